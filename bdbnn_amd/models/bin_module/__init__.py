from . import binarized_modules
