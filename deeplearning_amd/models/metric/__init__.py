from . import bdb  # noqa: F401
from .bdb import TripletLoss, hard_example_mining, pairwise_dist  # noqa: F401
from .evaluator import cmc_map  # noqa: F401
from .lovasz import lovasz_hinge, lovasz_softmax  # noqa: F401
