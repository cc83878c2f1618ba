from fengshen_amd.models.model_utils import (  # noqa: F401
    add_module_args,
    configure_optimizers,
    get_total_steps,
)
