from .rng import derive_seed, np_rng, torch_gen, sample_agents
from .logging import build_writer, print_exp_details, run_name
from .checkpoint import save_checkpoint, load_checkpoint
from .evaluation import get_loss_n_accuracy, materialize_eval_set

__all__ = ['derive_seed', 'np_rng', 'torch_gen', 'sample_agents',
           'build_writer', 'print_exp_details', 'run_name',
           'save_checkpoint', 'load_checkpoint', 'get_loss_n_accuracy',
           'materialize_eval_set']
