from .flash import flash_res_lse, hip_available, local_attention
from .reference import attention_reference

__all__ = ["flash_res_lse", "hip_available", "local_attention", "attention_reference"]
