from dlrover_amd.models.llama import LlamaConfig, LlamaForCausalLM  # noqa: F401
from dlrover_amd.models.nanogpt import GPTConfig, NanoGPT  # noqa: F401
