from .plots import ewma_vectorized, plot_run  # noqa: F401
