from .constants import *
