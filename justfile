# justfile — developer entry points (parity with the reference's justfile:
# run / querytest / test / test-all / docker builds; kind-e2e is replaced by
# the hermetic fake-apiserver e2e, which needs no cluster).

# dry-run the culler against a local/port-forwarded Prometheus, notebooks
# only, 48h idle window (mirrors the reference's `just run`)
run prometheus_url="http://localhost:9090":
    ./bin/gpu-pruner -e n -t 2880 --run-mode=dry-run --prometheus-url={{prometheus_url}}

# raw PromQL debug query
querytest query prometheus_url="http://localhost:9090":
    ./bin/querytest '{{query}}' {{prometheus_url}}

build:
    make -C native -j4 all

test: build
    make -C native unit
    python3 -m pytest tests -q -m "not gpu"

test-all: test
    make -C native asan
    make -C native tsan
    make -C native tsan-bin
    python3 -m pytest tests/test_tsan_binary.py -q

# GPU tier (requires an MI355X; the driver runs this via gpurun)
test-gpu:
    python3 -m pytest tests -q -m gpu

# real-apiserver tier: point KUBEBUILDER_ASSETS at a dir containing etcd +
# kube-apiserver (setup-envtest use -p path); skips cleanly otherwise
test-envtest:
    python3 -m pytest tests/test_envtest_e2e.py -q -rs

bench:
    python3 bench.py --steps 20 --warmup 3

build-docker:
    docker build -f deploy/Dockerfile -t gpu-pruner-amd:latest .
    docker build -f deploy/Dockerfile.exporter -t mi355-exporter:latest .

serve-exporter:
    ./bin/mi355-exporter --port 9400 --interval 1000
