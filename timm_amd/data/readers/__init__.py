from .reader_factory import create_reader
from .img_extensions import get_img_extensions, is_img_extension
