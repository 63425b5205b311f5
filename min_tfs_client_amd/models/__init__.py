"""Model families servable by the loopback PredictionService.

The reference serves TF SavedModels out of process; this framework serves
PyTorch-ROCm modules in process (random-init synthetic weights — no network
for checkpoints). Families match the BASELINE configs: ResNet-50 (config 2)
and BERT-base (config 3), plus the identity fixture model.
"""
from .resnet import resnet50, resnet50_servable  # noqa: F401
from .bert import bert_base, bert_servable  # noqa: F401
