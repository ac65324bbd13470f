# oracle/ — TEST INFRASTRUCTURE ONLY.
#
# CPU restatement of the reference hot-path arithmetic (Oobleck's fx-sharded
# GPT-2 layers: /root/reference/oobleck/module/sharding.py:12-47 layer
# boundaries, executed by /root/reference/oobleck/execution/pipeline.py:169-244).
#
# Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
# import this package, and only as the CHECKER / reported CPU baseline —
# never as the shipped compute path.  The product path (oobleck_amd) must
# fail loudly when the HIP extension is missing; it never falls back here.
from .gpt2_oracle import (  # noqa: F401
    OracleConfig,
    init_layer_params,
    layer_param_numel,
    layer_forward,
    model_forward,
    model_forward_backward,
    adamw_step,
)
