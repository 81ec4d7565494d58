"""Contract constants — parity with the reference's pkg/utils/consts
(SURVEY.md §8 ports/paths table)."""

PORT_INFERENCE = 5000            # OpenAI API (consts.go:140-141)
PORT_DECODE_ENGINE = 5001        # decode role engine behind router sidecar
PORT_KV_EVENTS = 5557            # KV event bus (consts.go:143-145)
PORT_RENDEZVOUS = 29500          # torchrun master (replaces Ray :6379)
PORT_TOKENIZER_SIDECAR = 8100
PORT_CONTROLLER_METRICS = 8080
PORT_CONTROLLER_PROBES = 8081

PATH_SHM = "/dev/shm"
PATH_CONFIG = "/mnt/config"
PATH_DATA = "/mnt/data"
PATH_ADAPTER = "/mnt/adapter"
PATH_WEIGHTS = "/workspace/weights"
NVME_STORAGE_CLASS = "kaito-local-nvme-disk"

ENV_INFERENCE_ROLE = "KAITO_INFERENCE_ROLE"
ENV_POD_INDEX = "POD_INDEX"
ENV_PROCESSOR = "KAITO_PROCESSOR"
ENV_ROCM_PATH = "ROCM_PATH"

BENCHMARK_CONFIG_TAG = "KAITO_BENCHMARK_CONFIG"
BENCHMARK_RESULT_TAG = "KAITO_BENCHMARK_RESULT"
