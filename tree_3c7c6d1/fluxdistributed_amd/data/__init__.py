from .loader import PrefetchLoader  # noqa: F401
from .synthetic import SyntheticBatcher  # noqa: F401
from .imagenet import minibatch, train_solutions, labels, makepaths  # noqa: F401
from .preprocess import preprocess, center_crop, resize_smallest_dimension  # noqa: F401
