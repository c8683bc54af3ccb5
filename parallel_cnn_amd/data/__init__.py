from .mnist import load_idx_images, load_idx_labels, load_mnist, synthetic_mnist  # noqa: F401
