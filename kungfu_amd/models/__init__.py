from kungfu_amd.models.slp import SLP  # noqa
from kungfu_amd.models.resnet import resnet50  # noqa
from kungfu_amd.models.bert import bert_base  # noqa
from kungfu_amd.models.vgg import vgg16  # noqa
from kungfu_amd.models.inception import inception_v3  # noqa
