from .bigfile_format import BigFile, FileMPI, Block, Dataset
