"""Pipeline runtime: a chained click group streaming task dicts through
generator operators.

Keeps the semantics of the reference's chunkflow/lib/flow.py: `main` is a
chain-mode click group (:44-59); each subcommand returns a function over the
task stream; the result callback threads one initial task through every
operator and drains the stream (:62-79); @operator/@generator mirror the
decorators at :82-105. A task is a plain dict carrying named chunks plus
'log': {'timer': {...}} and optionally 'bbox'/'bbox_index'.
"""
from functools import update_wrapper, wraps

import click

from .cartesian import Cartesian

# global pipeline state (reference lib/flow.py:20)
state = {'operators': {}}
DEFAULT_CHUNK_NAME = 'chunk'


def get_initial_task() -> dict:
    return {'log': {'timer': {}}}


def default_none(ctx, _, value):
    """click returns () for unset nargs=3 options; normalize to None
    (reference lib/flow.py:30-39)."""
    return value if value else None


class CartesianParamType(click.ParamType):
    name = 'Cartesian'

    def convert(self, value, param, ctx):
        assert len(value) == 3
        return Cartesian.from_collection(value)


CartesianParam = CartesianParamType()


@click.group(chain=True)
@click.option('--mip', '-m', type=click.INT, default=0,
              help='default mip level of chunks.')
@click.option('--dry-run/--real-run', default=False,
              help='dry run or real run. default is real run.')
@click.option('--verbose/--quiet', default=False,
              help='show more information or not. default is False.')
def main(mip, dry_run, verbose):
    """Compose operators and create your own pipeline."""
    state['mip'] = mip
    state['dry_run'] = dry_run
    state['verbose'] = verbose
    if dry_run:
        print('\nYou are using dry-run mode, will not do the work!')


@main.result_callback()
def process_commands(operators, mip, dry_run, verbose):
    stream = [get_initial_task()]
    for op in operators:
        stream = op(stream)
    for _ in stream:
        pass


def operator(func):
    """Subcommand body -> a function over the task stream."""
    @wraps(func)
    def wrapper(*args, **kwargs):
        def op(stream):
            return func(stream, *args, **kwargs)
        return op
    return wrapper


def generator(func):
    """Like operator, but the body yields fresh tasks and ignores the
    incoming stream contents."""
    @operator
    def new_func(stream, *args, **kwargs):
        for item in func(*args, **kwargs):
            yield item
    return update_wrapper(new_func, func)
