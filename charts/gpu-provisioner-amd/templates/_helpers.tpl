{{- define "gpu-provisioner-amd.name" -}}
{{- default .Chart.Name .Values.nameOverride | trunc 63 | trimSuffix "-" -}}
{{- end -}}

{{- define "gpu-provisioner-amd.fullname" -}}
{{- if .Values.fullnameOverride -}}
{{- .Values.fullnameOverride | trunc 63 | trimSuffix "-" -}}
{{- else -}}
{{- printf "%s" (include "gpu-provisioner-amd.name" .) -}}
{{- end -}}
{{- end -}}

{{- define "gpu-provisioner-amd.labels" -}}
helm.sh/chart: {{ printf "%s-%s" .Chart.Name .Chart.Version }}
app.kubernetes.io/name: {{ include "gpu-provisioner-amd.name" . }}
app.kubernetes.io/instance: {{ .Release.Name }}
app.kubernetes.io/version: {{ .Chart.AppVersion }}
app.kubernetes.io/managed-by: {{ .Release.Service }}
{{- end -}}

{{- define "gpu-provisioner-amd.selectorLabels" -}}
app.kubernetes.io/name: {{ include "gpu-provisioner-amd.name" . }}
app.kubernetes.io/instance: {{ .Release.Name }}
{{- end -}}

{{- define "gpu-provisioner-amd.serviceAccountName" -}}
{{- if .Values.serviceAccount.create -}}
{{- default (include "gpu-provisioner-amd.fullname" .) .Values.serviceAccount.name -}}
{{- else -}}
{{- default "default" .Values.serviceAccount.name -}}
{{- end -}}
{{- end -}}
