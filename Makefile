# gpu-provisioner-amd — target surface mirroring the reference Makefile
# (build, unit-test, e2etests, docker-build, az-* setup) adapted to the
# Python+HIP toolchain.

VERSION ?= 0.1.0
IMAGE ?= ghcr.io/kaito-project/gpu-provisioner-amd:$(VERSION)
PYTHON ?= python
HIPCC ?= hipcc
OFFLOAD_ARCH ?= gfx950

AZURE_SUBSCRIPTION_ID ?=
AZURE_RESOURCE_GROUP ?= gpu-provisioner-amd-rg
AZURE_CLUSTER_NAME ?= gpu-provisioner-amd-aks
AZURE_LOCATION ?= eastus2
IDENTITY_NAME ?= gpu-provisioner-amd-id

.PHONY: all build nodeagent unit-test gpu-test e2etests bench lint lint-full clean \
        docker-build docker-build-multiarch helm-template \
        az-mkrg az-mkaks az-identity az-federated-credential az-patch-helm

all: build

build: nodeagent

nodeagent: gpu_provisioner_amd/_native/libmi355x_nodeagent.so

gpu_provisioner_amd/_native/libmi355x_nodeagent.so: nodeagent/agent.hip
	mkdir -p gpu_provisioner_amd/_native
	$(HIPCC) --offload-arch=$(OFFLOAD_ARCH) -O3 -shared -fPIC $< -o $@

unit-test:
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

gpu-test: nodeagent
	$(PYTHON) -m pytest tests/ -q -m gpu

# the e2e suite: in-process backend, then the same 8 reference specs over
# the production HTTP transport (envtest-lite REST server)
e2etests:
	$(PYTHON) -m pytest tests/test_e2e_suite.py -q
	E2E_TRANSPORT=http $(PYTHON) -m pytest tests/test_e2e_suite.py -q

# live-cluster backend: requires E2E_LIVE=1 + KUBECONFIG (fails loudly without)
e2etests-live:
	E2E_LIVE=1 $(PYTHON) -m pytest tests/test_e2e_suite.py -q --timeout 3600

# reliability tiers: chaos (CHAOS_SEED=N sweeps interleavings), crash-restart,
# HA failover, workqueue/informer model checks
chaos-test:
	$(PYTHON) -m pytest tests/test_chaos.py tests/test_crash_restart.py \
	  tests/test_ha_failover.py tests/test_workqueue_model.py \
	  tests/test_informer_consistency.py -q

bench:
	$(PYTHON) bench.py --steps 10 --warmup 3

# offline tier: self-contained AST linter + bytecode compile; CI adds
# ruff + mypy on top (.github/workflows/lint.yml). `make lint-full` runs
# those too when they are installed locally.
lint:
	$(PYTHON) hack/lint.py
	$(PYTHON) -m compileall -q gpu_provisioner_amd tests hack bench.py

lint-full: lint
	@command -v ruff >/dev/null && ruff check . || echo "ruff not installed; skipped (CI runs it)"
	@command -v mypy >/dev/null && mypy gpu_provisioner_amd || echo "mypy not installed; skipped (CI runs it)"

docker-build:
	docker build --build-arg VERSION=$(VERSION) -t $(IMAGE) .

# multi-arch (reference Makefile:127-160 docker-buildx): PUSH=--push to publish
PLATFORMS ?= linux/amd64,linux/arm64
PUSH ?=
docker-build-multiarch:
	docker buildx build --platform $(PLATFORMS) $(PUSH) \
	  --build-arg VERSION=$(VERSION) -t $(IMAGE) .

helm-template:
	helm template gpu-provisioner-amd charts/gpu-provisioner-amd \
	  --namespace gpu-provisioner \
	  --set settings.azure.clusterName=$(AZURE_CLUSTER_NAME)

clean:
	rm -rf gpu_provisioner_amd/_native/*.so build/ .pytest_cache

# --- Azure cluster/identity setup (reference Makefile:62-122) ---------------

az-mkrg:
	az group create --name $(AZURE_RESOURCE_GROUP) --location $(AZURE_LOCATION)

az-mkaks:
	az aks create --name $(AZURE_CLUSTER_NAME) --resource-group $(AZURE_RESOURCE_GROUP) \
	  --node-count 1 --generate-ssh-keys --enable-oidc-issuer --enable-workload-identity

az-identity:
	az identity create --name $(IDENTITY_NAME) --resource-group $(AZURE_RESOURCE_GROUP)
	az role assignment create \
	  --assignee $$(az identity show --name $(IDENTITY_NAME) --resource-group $(AZURE_RESOURCE_GROUP) --query principalId -o tsv) \
	  --role "Contributor" \
	  --scope /subscriptions/$(AZURE_SUBSCRIPTION_ID)/resourceGroups/$(AZURE_RESOURCE_GROUP)

az-federated-credential:
	az identity federated-credential create --name gpu-provisioner-amd \
	  --identity-name $(IDENTITY_NAME) --resource-group $(AZURE_RESOURCE_GROUP) \
	  --issuer $$(az aks show --name $(AZURE_CLUSTER_NAME) --resource-group $(AZURE_RESOURCE_GROUP) --query oidcIssuerProfile.issuerUrl -o tsv) \
	  --subject system:serviceaccount:gpu-provisioner:gpu-provisioner-amd \
	  --audiences api://AzureADTokenExchange

az-patch-helm:
	./hack/deploy/configure-helm-values.sh $(AZURE_CLUSTER_NAME) $(AZURE_RESOURCE_GROUP) $(IDENTITY_NAME)
