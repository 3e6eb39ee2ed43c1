"""MAE self-supervised pretraining (ViT-B/16 encoder on visible patches,
8-block decoder, normalized-pixel MSE).  Reference capability: projects/MAE."""

from libai_amd.config import LazyCall
from libai_amd.data import build_image_train_loader
from libai_amd.data.datasets import SyntheticImageDataset
from libai_amd.models import MAEForPreTraining
from libai_amd.scheduler import WarmupCosineLR

from .common.optim import optim  # noqa: F401
from .common.train import train

model = LazyCall(MAEForPreTraining)(
    img_size=224,
    patch_size=16,
    embed_dim=768,
    depth=12,
    num_heads=12,
    decoder_embed_dim=512,
    decoder_depth=8,
    decoder_num_heads=16,
    mask_ratio=0.75,
    norm_pix_loss=True,
)

dataloader = dict(
    train=LazyCall(build_image_train_loader)(
        dataset=LazyCall(SyntheticImageDataset)(
            size=65536,
            img_size=224,
            num_classes=1000,
        ),
        train_batch_size=64,
        num_workers=4,
    ),
)

optim.lr = 1.5e-4
train.scheduler = LazyCall(WarmupCosineLR)(
    max_iter=2000,
    warmup_iter=200,
    warmup_factor=0.001,
    alpha=0.0,
)

train.update(
    output_dir="./output/mae_pretrain",
    train_micro_batch_size=64,
    train_iter=2000,
    log_period=10,
    amp=dict(enabled=True),
    evaluation=dict(enabled=False, eval_period=0),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
    ),
)
