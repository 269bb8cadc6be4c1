"""Plotting helpers for the examples (capability parity with reference
examples/plot.py: trimesh / vertex-scalar rendering and convergence-history
plots).  matplotlib only; safe to import headless (Agg backend).
"""
import math

import matplotlib

matplotlib.use("Agg")
import matplotlib.pyplot as plt  # noqa: E402
import numpy as np  # noqa: E402


def trimesh(vertices, indices, labels=False, ax=None):
    """Draw a 2-D triangle mesh; optionally label vertex ids."""
    from matplotlib import collections

    vertices = np.asarray(vertices)
    indices = np.asarray(indices)
    tris = vertices[indices.ravel(), :].reshape(
        (indices.shape[0], indices.shape[1], 2))
    col = collections.PolyCollection(
        tris, lw=1, edgecolor="black", facecolor="gray", alpha=0.5)
    if ax is None:
        _, ax = plt.subplots()
    ax.add_collection(col, autolim=True)
    ax.autoscale_view()
    if labels:
        for i, (x, y) in enumerate(vertices):
            ax.annotate(str(i), (x, y))
    return ax


def vertex_scalar(vertices, values, ax=None, cmap="viridis", s=4):
    """Scatter a per-vertex scalar field over 2-D vertex positions."""
    vertices = np.asarray(vertices)
    values = np.asarray(values)
    if ax is None:
        _, ax = plt.subplots()
    sc = ax.scatter(vertices[:, 0], vertices[:, 1], c=values, cmap=cmap, s=s)
    plt.colorbar(sc, ax=ax)
    return ax


def grid_scalar(values, nx=None, ax=None, cmap="viridis"):
    """Render a flattened nx*ny grid field (e.g. a Poisson solution)."""
    values = np.asarray(values)
    if nx is None:
        nx = int(round(math.sqrt(values.size)))
    img = values.reshape(-1, nx)
    if ax is None:
        _, ax = plt.subplots()
    im = ax.imshow(img, origin="lower", cmap=cmap)
    plt.colorbar(im, ax=ax)
    return ax


def convergence_history(residuals, labels=None, ax=None, fname=None):
    """Semilog plot of one or more residual histories."""
    if ax is None:
        _, ax = plt.subplots()
    rs = residuals if isinstance(residuals, (list, tuple)) and \
        hasattr(residuals[0], "__len__") else [residuals]
    for i, r in enumerate(rs):
        lbl = labels[i] if labels else f"run {i}"
        ax.semilogy(np.asarray(r), label=lbl)
    ax.set_xlabel("iteration")
    ax.set_ylabel("residual norm")
    ax.legend()
    if fname:
        ax.figure.savefig(fname, bbox_inches="tight", dpi=120)
    return ax
