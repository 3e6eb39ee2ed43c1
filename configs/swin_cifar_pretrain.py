"""Swin-T image classification on synthetic images.

Reference recipe: configs/swin_cifar100.py (SwinTransformer, mixup, AdamW).
"""

from libai_amd.config import LazyCall
from libai_amd.data import build_image_train_loader
from libai_amd.data.datasets import SyntheticImageDataset
from libai_amd.data.mixup import Mixup
from libai_amd.models import SwinTransformer

from .common.optim import optim
from .common.train import train

model = LazyCall(SwinTransformer)(
    img_size=224,
    patch_size=4,
    embed_dim=96,
    depths=(2, 2, 6, 2),
    num_heads=(3, 6, 12, 24),
    window_size=7,
    num_classes=100,
    drop_path_rate=0.2,
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
    output_dir="./output/swin_pretrain",
    train_micro_batch_size=64,
    train_iter=1000,
    log_period=10,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=12,
    ),
)
