"""Real-data acceptance: meta-train MAML++ on actual Omniglot episodes
(compact 28x28 npz shipped in-repo) on the GPU and check few-shot val
accuracy.  This is the accuracy-parity smoke for the full stack: real
episode sampler + bf16 HIP kernels + second-order + MSL + LSLR."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from howtotrainyourmamlpytorch_amd.config import get_args
from howtotrainyourmamlpytorch_amd.data import MetaLearningSystemDataLoader
from howtotrainyourmamlpytorch_amd.meta.engine import MAMLFewShotClassifier


def test_omniglot_5way_1shot_quick_training():
    args = get_args([
        "--dataset_name", "omniglot_dataset",
        "--dataset_path", "datasets/omniglot_28x28.npz",
        "--batch_size", "16",
        "--num_classes_per_set", "5",
        "--num_samples_per_class", "1",
        "--num_target_samples", "1",
        "--image_height", "28", "--image_width", "28", "--image_channels", "1",
        "--cnn_num_filters", "64",
        "--number_of_training_steps_per_iter", "5",
        "--number_of_evaluation_steps_per_iter", "5",
        "--multi_step_loss_num_epochs", "10",
        "--total_epochs", "100", "--total_iter_per_epoch", "500",
        "--meta_learning_rate", "0.001",
        "--num_dataprovider_workers", "2",
        "--seed", "104",
    ])
    device = torch.device("cuda", 0)
    model = MAMLFewShotClassifier(im_shape=(2, 1, 28, 28), device=device, args=args)
    loader = MetaLearningSystemDataLoader(args)

    accs = []
    for i, batch in enumerate(loader.get_train_batches(300, augment_images=True)):
        losses, _ = model.run_train_iter(batch, epoch=0)
        accs.append(losses["accuracy"])
    early = sum(accs[:20]) / 20
    late = sum(accs[-20:]) / 20

    val_accs = []
    for batch in loader.get_val_batches(8):
        vl, _ = model.run_validation_iter(batch)
        val_accs.append(vl["accuracy"])
    val_acc = sum(val_accs) / len(val_accs)
    print(f"omniglot 5w1s: train early={early:.3f} late={late:.3f} val={val_acc:.3f}")
    # 5-way chance = 0.2; 300 iters of MAML++ on Omniglot must be well
    # into learning (full runs reach ~99%)
    assert late > 0.75, f"train acc too low: early={early:.3f} late={late:.3f}"
    assert val_acc > 0.70, f"val acc too low: {val_acc:.3f}"
