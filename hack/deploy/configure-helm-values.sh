#!/usr/bin/env bash
# Render gpu-provisioner-amd-values.yaml from the live AKS cluster + managed
# identity (reference hack/deploy/configure-helm-values.sh:1-35).
set -euo pipefail

CLUSTER_NAME=${1:?usage: $0 <cluster-name> <resource-group> <identity-name>}
RESOURCE_GROUP=${2:?usage: $0 <cluster-name> <resource-group> <identity-name>}
IDENTITY_NAME=${3:?usage: $0 <cluster-name> <resource-group> <identity-name>}

SUBSCRIPTION_ID=$(az account show --query id -o tsv)
TENANT_ID=$(az account show --query tenantId -o tsv)
LOCATION=$(az aks show --name "$CLUSTER_NAME" --resource-group "$RESOURCE_GROUP" --query location -o tsv)
IDENTITY_CLIENT_ID=$(az identity show --name "$IDENTITY_NAME" --resource-group "$RESOURCE_GROUP" --query clientId -o tsv)

cat > gpu-provisioner-amd-values.yaml <<EOF
settings:
  azure:
    location: "${LOCATION}"
    resourceGroup: "${RESOURCE_GROUP}"
    subscriptionID: "${SUBSCRIPTION_ID}"
    tenantID: "${TENANT_ID}"
    clusterName: "${CLUSTER_NAME}"
    clientID: "${IDENTITY_CLIENT_ID}"
  deploymentMode: self-hosted
serviceAccount:
  annotations:
    azure.workload.identity/client-id: "${IDENTITY_CLIENT_ID}"
    azure.workload.identity/tenant-id: "${TENANT_ID}"
EOF

echo "wrote gpu-provisioner-amd-values.yaml"
