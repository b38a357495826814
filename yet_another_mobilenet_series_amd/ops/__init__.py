from .backend import (  # noqa: F401
    engine_mode,
    extension_available,
    load_extension,
    set_engine_mode,
    use_hip,
)
from .functional import (  # noqa: F401
    ACT_HSWISH,
    ACT_IDENTITY,
    ACT_RELU,
    ACT_RELU6,
    conv_bn_act,
    global_avg_pool,
    label_smooth_ce,
    linear,
    squeeze_excite,
)
