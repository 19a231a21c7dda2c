"""Drop-in import shim for the reference's utils package."""
from . import utils
from . import KD_loss
