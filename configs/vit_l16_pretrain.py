"""ViT-L/16 ImageNet-shape synthetic (BASELINE config #5: TP=2 PP=4)."""

from libai_amd.config import ConfigDict, LazyCall
from libai_amd.data import build_image_train_loader
from libai_amd.data.datasets import SyntheticImageDataset
from libai_amd.models import VisionTransformer

from .common.optim import optim
from .common.train import train

vit_cfg = ConfigDict(
    img_size=224,
    patch_size=16,
    embed_dim=1024,
    depth=24,
    num_heads=16,
    mlp_ratio=4.0,
    num_classes=1000,
)

model = LazyCall(VisionTransformer)(cfg=vit_cfg)

dataloader = dict(
    train=LazyCall(build_image_train_loader)(
        dataset=LazyCall(SyntheticImageDataset)(img_size=224, num_classes=1000,
                                                size=65536),
        train_batch_size=32,
        num_workers=2,
    ),
)

train.update(
    output_dir="./output/vit_l16_pretrain",
    train_micro_batch_size=32,
    num_accumulation_steps=4,
    train_iter=1000,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=1,
        tensor_parallel_size=2,
        pipeline_parallel_size=4,
        pipeline_num_layers=vit_cfg.depth,
    ),
)
