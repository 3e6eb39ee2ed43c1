"""ConvNeXt-T image classification on synthetic images.

Reference recipe: projects/ConvNeXT (ConvNeXt on the library; DP training,
mixup, AdamW — same training harness as the Swin recipe).
"""

from libai_amd.config import LazyCall
from libai_amd.data import build_image_train_loader
from libai_amd.data.datasets import SyntheticImageDataset
from libai_amd.data.mixup import Mixup
from libai_amd.models import ConvNeXt

from .common.optim import optim
from .common.train import train

model = LazyCall(ConvNeXt)(
    img_size=224,
    num_classes=100,
    depths=(3, 3, 9, 3),
    dims=(96, 192, 384, 768),
    drop_path_rate=0.1,
)

dataloader = dict(
    train=LazyCall(build_image_train_loader)(
        dataset=LazyCall(SyntheticImageDataset)(size=8192, img_size=224,
                                                num_classes=100),
        train_batch_size=64,
        num_workers=2,
        mixup_func=LazyCall(Mixup)(mixup_alpha=0.8, cutmix_alpha=1.0,
                                   label_smoothing=0.1, num_classes=100),
    ),
)

train.update(
    output_dir="./output/convnext_pretrain",
    train_micro_batch_size=64,
    train_iter=1000,
    log_period=10,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=18,
    ),
)
