from .fftpower import FFTPower, FFTBase, project_to_basis
