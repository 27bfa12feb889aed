from .gpt2 import GPT2Config, GPT2ForCausalLM
from .llama import LlamaConfig, LlamaForCausalLM
from .lora import (FrozenLinear, LoRALinearModule, load_adapter,
                   lora_state_dict, save_adapter)

MODEL_REGISTRY = {
    "llama2-7b": lambda **kw: LlamaForCausalLM(LlamaConfig.llama2_7b(), **kw),
    "llama2-13b": lambda **kw: LlamaForCausalLM(LlamaConfig.llama2_13b(), **kw),
    "gpt2-small": lambda **kw: GPT2ForCausalLM(GPT2Config.small(), **kw),
}
