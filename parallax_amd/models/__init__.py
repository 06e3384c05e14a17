from parallax_amd.models.simple import build_simple_graph
from parallax_amd.models.lm1b import LM1BModel, build_lm1b_graph
from parallax_amd.models.resnet import ResNet, build_resnet_graph
from parallax_amd.models.cnn import MODEL_REGISTRY, build_cnn_graph
from parallax_amd.models.nmt import (NMTTransformer, GNMTAttention,
                                     build_nmt_graph)
from parallax_amd.models.skip_thoughts import (SkipThoughts,
                                               build_skip_thoughts_graph)
