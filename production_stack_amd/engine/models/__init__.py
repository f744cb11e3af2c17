from production_stack_amd.engine.models.llama import LlamaForCausalLM

__all__ = ["LlamaForCausalLM"]
