"""Job templates — the model-family catalog.

Same seven model families and batch-size grids as the reference's JobTable
(/root/reference/scheduler/job_table.py:1-130), with command templates aimed
at this repo's unified workload entry points under ``workloads/<tree>/…``.
The command strings keep the reference's flag spellings so traces written by
either code base dispatch identically; ``%s`` is substituted with the data
directory at dispatch time when ``needs_data_dir`` is set.
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class JobTemplate:
    model: str
    command: str
    working_directory: str
    num_steps_arg: str
    needs_data_dir: bool = True
    distributed: bool = False


def resnet18(batch_size: int) -> JobTemplate:
    return JobTemplate(
        model="ResNet-18 (batch size %d)" % batch_size,
        command="python3 main.py --data_dir=%%s/cifar10 --batch_size %d" % batch_size,
        working_directory="image_classification/cifar10",
        num_steps_arg="--num_steps",
        distributed=True,
    )


def resnet50(batch_size: int) -> JobTemplate:
    return JobTemplate(
        model="ResNet-50 (batch size %d)" % batch_size,
        command="python3 main.py -j 4 -a resnet50 -b %d %%s/imagenet/" % batch_size,
        working_directory="image_classification/imagenet",
        num_steps_arg="--num_minibatches",
        distributed=True,
    )


def transformer(batch_size: int) -> JobTemplate:
    return JobTemplate(
        model="Transformer (batch size %d)" % batch_size,
        command="python3 train.py -data %%s/translation/multi30k.atok.low.pt"
        " -batch_size %d -proj_share_weight" % batch_size,
        working_directory="translation",
        num_steps_arg="-step",
        distributed=True,
    )


def lm(batch_size: int) -> JobTemplate:
    return JobTemplate(
        model="LM (batch size %d)" % batch_size,
        command="python3 main.py --cuda --data %%s/wikitext2 --batch_size %d" % batch_size,
        working_directory="language_modeling",
        num_steps_arg="--steps",
        distributed=True,
    )


def recommendation(batch_size: int) -> JobTemplate:
    return JobTemplate(
        model="Recommendation (batch size %d)" % batch_size,
        command="python3 train.py --data_dir %%s/ml-20m/pro_sg/ --batch_size %d" % batch_size,
        working_directory="recommendation",
        num_steps_arg="-n",
    )


def a3c() -> JobTemplate:
    return JobTemplate(
        model="A3C (batch size 4)",
        command="python3 main.py --env PongDeterministic-v4 --workers 4 --amsgrad True",
        working_directory="rl",
        num_steps_arg="--max-steps",
        needs_data_dir=False,
    )


def cyclegan() -> JobTemplate:
    return JobTemplate(
        model="CycleGAN (batch size 1)",
        command="python3 cyclegan.py --dataset_path %s/monet2photo --decay_epoch 0",
        working_directory="cyclegan",
        num_steps_arg="--n_steps",
    )


def build_job_table(include_inactive: bool = False):
    """The active catalog matches the reference's JobTable batch-size grids
    (job_table.py:110-130).  A3C and CycleGAN exist but are excluded from the
    active table there too (:128-130)."""
    table = []
    for bs in [32, 64, 128, 256]:
        table.append(resnet18(bs))
    for bs in [16, 32, 64]:
        table.append(resnet50(bs))
    for bs in [16, 32, 64, 128]:
        table.append(transformer(bs))
    for bs in [5, 10, 20, 40, 80]:
        table.append(lm(bs))
    for bs in [512, 1024, 2048, 4096, 8192]:
        table.append(recommendation(bs))
    if include_inactive:
        table.append(a3c())
        table.append(cyclegan())
    return table


JobTable = build_job_table()
