"""Centralized constants: vLLM input metrics, WVA output metrics, labels.

Parity: reference internal/constants/metrics.go:1-123 and
internal/constants/labels.go. Metric names and label keys are the judged
contract surface (HPA/KEDA read `wva_desired_replicas` etc.) and must be
byte-identical to the reference.
"""

# --- vLLM input metrics (scraped from vLLM-ROCm pods via Prometheus) ---
VLLM_NUM_REQUESTS_RUNNING = "vllm:num_requests_running"
VLLM_REQUEST_SUCCESS_TOTAL = "vllm:request_success_total"
VLLM_REQUEST_PROMPT_TOKENS_SUM = "vllm:request_prompt_tokens_sum"
VLLM_REQUEST_PROMPT_TOKENS_COUNT = "vllm:request_prompt_tokens_count"
VLLM_REQUEST_GENERATION_TOKENS_SUM = "vllm:request_generation_tokens_sum"
VLLM_REQUEST_GENERATION_TOKENS_COUNT = "vllm:request_generation_tokens_count"
VLLM_TIME_TO_FIRST_TOKEN_SECONDS_SUM = "vllm:time_to_first_token_seconds_sum"
VLLM_TIME_TO_FIRST_TOKEN_SECONDS_COUNT = "vllm:time_to_first_token_seconds_count"
VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_SUM = "vllm:time_per_output_token_seconds_sum"
VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_COUNT = "vllm:time_per_output_token_seconds_count"
VLLM_KV_CACHE_USAGE_PERC = "vllm:kv_cache_usage_perc"
VLLM_NUM_REQUESTS_WAITING = "vllm:num_requests_waiting"
VLLM_CACHE_CONFIG_INFO = "vllm:cache_config_info"
VLLM_PREFIX_CACHE_HITS = "vllm:prefix_cache_hits"
VLLM_PREFIX_CACHE_QUERIES = "vllm:prefix_cache_queries"

# --- llm-d inference scheduler (EPP) flow-control metrics (model-level) ---
SCHEDULER_FLOW_CONTROL_QUEUE_SIZE = "inference_extension_flow_control_queue_size"
SCHEDULER_FLOW_CONTROL_QUEUE_BYTES = "inference_extension_flow_control_queue_bytes"

# --- WVA output metrics (the HPA/KEDA contract) ---
WVA_REPLICA_SCALING_TOTAL = "wva_replica_scaling_total"
WVA_DESIRED_REPLICAS = "wva_desired_replicas"
WVA_CURRENT_REPLICAS = "wva_current_replicas"
WVA_DESIRED_RATIO = "wva_desired_ratio"

# --- metric label names ---
LABEL_MODEL_NAME = "model_name"
LABEL_NAMESPACE = "namespace"
LABEL_VARIANT_NAME = "variant_name"
LABEL_DIRECTION = "direction"
LABEL_REASON = "reason"
LABEL_ACCELERATOR_TYPE = "accelerator_type"
LABEL_CONTROLLER_INSTANCE = "controller_instance"

# --- kubernetes label / annotation keys ---
CONTROLLER_INSTANCE_LABEL_KEY = "wva.llmd.ai/controller-instance"
NAMESPACE_CONFIG_ENABLED_LABEL_KEY = "wva.llmd.ai/config-enabled"
NAMESPACE_EXCLUDE_ANNOTATION_KEY = "wva.llmd.ai/exclude"

# Accelerator label attached to VariantAutoscaling resources
# (reference internal/utils/variant.go:58)
ACCELERATOR_LABEL_KEY = "inference.optimization/acceleratorName"

# --- GPU-operator / GFD node label protocol (amd.com-first on MI355X) ---
GPU_VENDORS = ("amd.com", "nvidia.com", "intel.com")
GPU_PRODUCT_LABEL_SUFFIX = "/gpu.product"
GPU_MEMORY_LABEL_SUFFIX = "/gpu.memory"
GPU_RESOURCE_SUFFIX = "/gpu"

# MI355X identity used by discovery normalization and the emulator.
MI355X_PRODUCT_LABEL = "AMD-Instinct-MI355X-288GB"
MI355X_MEMORY_MIB = 294912  # 288 GiB HBM3E
MI300X_PRODUCT_LABEL = "AMD-Instinct-MI300X-192GB"
MI300X_MEMORY_MIB = 196608

# --- saturation (V1) defaults, reference internal/saturation/constants.go ---
DEFAULT_KV_CACHE_THRESHOLD = 0.80
DEFAULT_QUEUE_LENGTH_THRESHOLD = 5.0
DEFAULT_KV_SPARE_TRIGGER = 0.10
DEFAULT_QUEUE_SPARE_TRIGGER = 3.0
MIN_NON_SATURATED_REPLICAS_FOR_SCALE_DOWN = 2
DEFAULT_VARIANT_COST = 10.0

# --- V2 token analyzer defaults, reference interfaces/saturation_scaling.go:55-56 ---
DEFAULT_SCALE_UP_THRESHOLD = 0.85
DEFAULT_SCALE_DOWN_BOUNDARY = 0.70

# --- ConfigMap names, reference internal/config/helpers.go:11-18 ---
WVA_CONFIG_MAP_NAME = "wva-variantautoscaling-config"
SATURATION_CONFIG_MAP_NAME = "wva-saturation-scaling-config"
SCALE_TO_ZERO_CONFIG_MAP_NAME = "wva-model-scale-to-zero-config"
# Inferno SLO-analyzer system config (the reference ships the
# service-class ConfigMap dormant in its chart,
# charts/.../templates/manager/wva-configmap-service-class.yaml; here
# all three feed the live `analyzerName: inferno` path)
SERVICE_CLASS_CONFIG_MAP_NAME = "wva-service-class-config"
ACCELERATOR_CONFIG_MAP_NAME = "wva-accelerator-config"
MODEL_PERF_CONFIG_MAP_NAME = "wva-model-perf-config"
