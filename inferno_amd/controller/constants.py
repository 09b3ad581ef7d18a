"""Metric-name constants — byte-identical to the reference
(internal/constants/metrics.go:7-79): the vLLM input series queried from
Prometheus and the inferno_* output series consumed by HPA/KEDA."""

# vLLM input metrics
VLLM_NUM_REQUEST_RUNNING = "vllm:num_requests_running"
VLLM_REQUEST_SUCCESS_TOTAL = "vllm:request_success_total"
VLLM_REQUEST_PROMPT_TOKENS_SUM = "vllm:request_prompt_tokens_sum"
VLLM_REQUEST_PROMPT_TOKENS_COUNT = "vllm:request_prompt_tokens_count"
VLLM_REQUEST_GENERATION_TOKENS_SUM = "vllm:request_generation_tokens_sum"
VLLM_REQUEST_GENERATION_TOKENS_COUNT = "vllm:request_generation_tokens_count"
VLLM_TIME_TO_FIRST_TOKEN_SECONDS_SUM = "vllm:time_to_first_token_seconds_sum"
VLLM_TIME_TO_FIRST_TOKEN_SECONDS_COUNT = "vllm:time_to_first_token_seconds_count"
VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_SUM = "vllm:time_per_output_token_seconds_sum"
VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_COUNT = "vllm:time_per_output_token_seconds_count"

# Inferno output metrics
INFERNO_REPLICA_SCALING_TOTAL = "inferno_replica_scaling_total"
INFERNO_DESIRED_REPLICAS = "inferno_desired_replicas"
INFERNO_CURRENT_REPLICAS = "inferno_current_replicas"
INFERNO_DESIRED_RATIO = "inferno_desired_ratio"

# label names
LABEL_MODEL_NAME = "model_name"
LABEL_NAMESPACE = "namespace"
LABEL_VARIANT_NAME = "variant_name"
LABEL_DIRECTION = "direction"
LABEL_REASON = "reason"
LABEL_ACCELERATOR_TYPE = "accelerator_type"
