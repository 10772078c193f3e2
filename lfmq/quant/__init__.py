from .rtn import RTNQuantizer
from .gptq import GPTQQuantizer, gptq_quantize_layer
from .awq import AWQQuantizer, search_scale, apply_scale
from .calib import CalibStream, InputCatcher
