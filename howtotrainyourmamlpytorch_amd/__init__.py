"""MI355X-native MAML / MAML++ few-shot meta-learning framework.

A from-scratch re-design of the capabilities of
AntreasAntoniou/HowToTrainYourMAMLPytorch (see /root/reference and SURVEY.md)
for AMD Instinct MI355X (CDNA4 / gfx950):

* the whole meta-batch of inner loops runs **task-batched** in one autograd
  graph over a flat fast-weight arena ``[tasks, P]`` (the reference loops
  tasks serially in Python, ``few_shot_learning_system.py:193``);
* the hot ops are hand-written HIP/CDNA4 kernels (MFMA 3x3 conv, fused
  per-step BN + leaky-ReLU, fused LSLR arena update, fused softmax-CE,
  fused Adam) with a pure-PyTorch reference path used on CPU and as the
  numerics oracle;
* multi-GPU runs are one process per GPU with a single flat RCCL
  all-reduce of meta-gradients over xGMI (the reference uses
  ``nn.DataParallel`` over the within-task image batch,
  ``few_shot_learning_system.py:77``).

The public CLI (``train_maml_system.py``), the ``experiment_config`` JSON
schema and the checkpoint layout stay compatible with the reference.
"""

__version__ = "0.1.0"
