from parallax_amd.models.simple import build_simple_graph
from parallax_amd.models.lm1b import LM1BModel, build_lm1b_graph
from parallax_amd.models.resnet import ResNet, build_resnet_graph
