from .qlinear import QLinear, Int8PerTensorLinear, ColumnParallelQLinear, RowParallelQLinear
from .llama import LlamaConfig, LlamaForCausalLM
from .opt import OPTConfig, OPTForCausalLM
from .kvcache import KVCache
