"""Main entrypoint for reproduction runs — phase CLI compatible with the
reference's reproduction.py (same --phase values, same case-study names,
same run-id semantics [-1 = all, 0..99]), plus non-interactive flags so it
can run headless (--case-study / --run / --yes / --eval-type).

Reference: /root/reference/reproduction.py:12-204.
"""

import logging
import os
from enum import Enum
from typing import Optional

import click
import typer

from simple_tip_amd.config import MAX_NUM_MODELS, OUTPUT_FOLDER


class ReproductionType(str, Enum):
    TRAINING = "training"
    TEST_PRIO = "test_prio"
    ACTIVE_LEARNING = "active_learning"
    EVAL = "evaluation"
    ACTIVATION_COLLECTION = "at_collection"


class CaseStudyType(str, Enum):
    MNIST = "mnist"
    CIFAR10 = "cifar10"
    FASHION_MNIST = "fmnist"
    IMDB = "imdb"
    CIFAR10_RESNET = "cifar10_resnet"


class EvalType(str, Enum):
    TEST_PRIO = "test_prio"
    ACTIVE_LEARNING = "active_learning"
    APFD_STATS = "test_prio_statistics"
    ACTIVE_STATS = "active_learning_statistics"


app = typer.Typer(add_completion=False)


def _run_eval(eval_type: str):
    from simple_tip_amd.results import apfd_table, active_table, correlation

    if eval_type == EvalType.TEST_PRIO.value:
        apfd_table.run()
    elif eval_type == EvalType.ACTIVE_LEARNING.value:
        active_table.run()
    elif eval_type == EvalType.APFD_STATS.value:
        correlation.run_apfd()
    elif eval_type == EvalType.ACTIVE_STATS.value:
        correlation.run_active()
    else:
        raise ValueError(f"Unknown eval type: {eval_type}")
    typer.echo(f"Done. Results under {OUTPUT_FOLDER}/results/")


def main(
    phase: ReproductionType = typer.Option(
        "evaluation", prompt="Please select the type of work to reproduce"
    ),
    case_study: Optional[CaseStudyType] = typer.Option(None),
    run: Optional[int] = typer.Option(
        None, help="Run id to reproduce; -1 for all runs [0..99]"
    ),
    eval_type: Optional[EvalType] = typer.Option(None),
    num_processes: int = typer.Option(0, help="0 = run inline"),
    yes: bool = typer.Option(False, "--yes", "-y", help="skip confirmations"),
):
    """Reproduce training / test-prio / active-learning / evaluation phases."""
    logging.basicConfig(level=logging.INFO)
    os.makedirs(OUTPUT_FOLDER, exist_ok=True)

    if phase == ReproductionType.EVAL:
        if eval_type is None:
            eval_type = typer.prompt(
                "Which outcome do you want to reproduce?",
                type=click.Choice([c.value for c in EvalType], case_sensitive=False),
            )
        _run_eval(eval_type.value if isinstance(eval_type, EvalType) else eval_type)
        return

    if case_study is None:
        case_study = typer.prompt(
            "Please enter the case study you want to run",
            type=click.Choice([c.value for c in CaseStudyType], case_sensitive=False),
        )
    cs_name = case_study.value if isinstance(case_study, CaseStudyType) else case_study
    if run is None:
        run = typer.prompt(
            "Please enter the run(s) you want to reproduce (-1 for all) [-1, 0-99]",
            type=int,
        )
    if run == -1:
        if not yes:
            typer.confirm(
                f"Reproduce all {MAX_NUM_MODELS} runs for {cs_name}? This may "
                "take a long time.",
                default=False,
                abort=True,
            )
        run_ids = list(range(MAX_NUM_MODELS))
    else:
        run_ids = [run]

    from simple_tip_amd.studies import get_case_study

    cs_runner = get_case_study(cs_name)
    if phase == ReproductionType.TRAINING:
        cs_runner.train(run_ids, num_processes=num_processes)
    elif phase == ReproductionType.TEST_PRIO:
        cs_runner.run_prio_eval(run_ids, num_processes=num_processes)
    elif phase == ReproductionType.ACTIVE_LEARNING:
        cs_runner.run_active_learning_eval(run_ids, num_processes=num_processes)
    elif phase == ReproductionType.ACTIVATION_COLLECTION:
        cs_runner.collect_activations(run_ids, num_processes=num_processes)
    else:
        raise ValueError(f"Unknown phase: {phase}")
    typer.echo("Done.")


if __name__ == "__main__":
    typer.run(main)
