from horovod_amd.models.resnet import (ResNet, resnet50, resnet101,  # noqa: F401
                                       resnet152)
from horovod_amd.models.mlp import MNISTNet  # noqa: F401
from horovod_amd.models.bert import (BertForPretraining, bert_base,  # noqa: F401
                                     bert_large)
