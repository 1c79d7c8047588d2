from .builders import (build_half_plus_two, build_mlp, build_resnet50,
                       build_bert, write_model_repo)  # noqa: F401
