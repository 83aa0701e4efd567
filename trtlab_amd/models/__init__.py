"""trtlab_amd.models — model-family IR builders (random-init weights,
synthetic-data protocol, matching the reference's models/README.md:4-8
'random weights ... only good for synthetic tests')."""
from trtlab_amd.models.bert import build_bert  # noqa: F401
from trtlab_amd.models.gpt2 import build_gpt2  # noqa: F401
from trtlab_amd.models.resnet import build_resnet  # noqa: F401
from trtlab_amd.models.llama import build_llama  # noqa: F401
from trtlab_amd.models.vit import build_vit  # noqa: F401
from trtlab_amd.models.densenet import build_densenet  # noqa: F401
from trtlab_amd.models.checkpoint import (  # noqa: F401
    build_gpt2_from_safetensors, build_llama_from_safetensors)
