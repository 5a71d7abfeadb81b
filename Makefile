# Dev workflow — the reference's Makefile roles (/root/reference/Makefile,
# /root/reference/acp/Makefile) for the MI355X-native build.

PY ?= python3
KIND_CLUSTER ?= acp-amd

.PHONY: help build test test-gpu bench serve fmt crds release-manifests \
        docker-build kind-up kind-down deploy-local-kind observability-up

help: ## list targets
	@grep -E '^[a-zA-Z_-]+:.*## ' $(MAKEFILE_LIST) | awk -F':.*## ' '{printf "  %-22s %s\n", $$1, $$2}'

build: ## compile the gfx950 HIP extension in-tree (cross-compiles without a GPU)
	$(PY) build_ext.py

test: ## CPU test suite (the driver's round check)
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu: ## kernel/engine tests vs fp32 oracles (needs an MI355X)
	$(PY) -m pytest tests -q -m gpu

bench: ## flagship agent-loop benchmark (BASELINE.json config 3)
	$(PY) bench.py --steps 10 --warmup 2

serve: ## control plane + engine + REST on :8082
	$(PY) -m agentcontrolplane_amd serve --auto-approve

crds: ## regenerate config/crd/bases + the release CRD manifest
	$(PY) tools/gen_crds.py

release-manifests: crds ## rebuild config/release/latest.yaml from parts
	$(PY) -c "parts=[open(p).read().strip() for p in ['config/release/latest-crd.yaml','config/rbac/service_account.yaml','config/rbac/role.yaml','config/rbac/role_binding.yaml','config/rbac/leader_election_role.yaml','config/rbac/leader_election_role_binding.yaml','config/manager/manager.yaml']]; open('config/release/latest.yaml','w').write('\n'.join(parts)+'\n')"

docker-build: ## controller image (ROCm base; bundles python for stdio MCP)
	docker build -t acp-amd-controller:latest -f deploy/Dockerfile .

kind-up: ## create a kind cluster (reference Makefile:44-120 flow)
	kind create cluster --name $(KIND_CLUSTER)

kind-down:
	kind delete cluster --name $(KIND_CLUSTER)

deploy-local-kind: docker-build ## build + load + install into kind
	kind load docker-image acp-amd-controller:latest --name $(KIND_CLUSTER)
	kubectl apply -k config/default

observability-up: ## collector -> Tempo + Prometheus + Grafana (acp-example role)
	docker compose -f observability/docker-compose.yaml up -d
