"""Visualization suite.

Replicates reference mrcnn/visualize.py (P18, ~500 LoC): display_images,
random_colors, apply_mask, display_instances, display_differences,
draw_rois, draw_box, display_top_masks, plot_precision_recall,
plot_overlaps, draw_boxes, display_table, display_weight_stats.
matplotlib-based; contour drawing uses a marching-squares implementation
in-repo (skimage is unavailable).
"""

from __future__ import annotations

import colorsys
import random

import numpy as np

from .utils.metrics import compute_matches


def _plt():
    import matplotlib
    matplotlib.use("Agg", force=False)
    import matplotlib.pyplot as plt
    return plt


def find_contours(mask: np.ndarray, level: float = 0.5):
    """Minimal marching-squares contour extraction (skimage stand-in).

    Returns a list of [N, (y, x)] arrays tracing boundaries of mask>level.
    Good enough for plotting; not topology-exact for all corner cases.
    """
    m = (mask > level).astype(np.uint8)
    H, W = m.shape
    segments = []
    # Boundary edges between binary cells.
    for y in range(H):
        for x in range(W):
            if not m[y, x]:
                continue
            if y == 0 or not m[y - 1, x]:
                segments.append(((y - 0.5, x - 0.5), (y - 0.5, x + 0.5)))
            if y == H - 1 or not m[y + 1, x]:
                segments.append(((y + 0.5, x - 0.5), (y + 0.5, x + 0.5)))
            if x == 0 or not m[y, x - 1]:
                segments.append(((y - 0.5, x - 0.5), (y + 0.5, x - 0.5)))
            if x == W - 1 or not m[y, x + 1]:
                segments.append(((y - 0.5, x + 0.5), (y + 0.5, x + 0.5)))
    if not segments:
        return []
    # Chain segments into polylines.
    from collections import defaultdict
    adj = defaultdict(list)
    for a, b in segments:
        adj[a].append(b)
        adj[b].append(a)
    visited = set()
    contours = []
    for a, b in segments:
        if (a, b) in visited or (b, a) in visited:
            continue
        path = [a, b]
        visited.add((a, b))
        cur, prev = b, a
        while True:
            nxts = [p for p in adj[cur]
                    if p != prev and (cur, p) not in visited and
                    (p, cur) not in visited]
            if not nxts:
                break
            nxt = nxts[0]
            visited.add((cur, nxt))
            path.append(nxt)
            prev, cur = cur, nxt
            if nxt == path[0]:
                break
        contours.append(np.array(path, dtype=np.float32))
    return contours


def random_colors(N, bright=True):
    """HSV-spaced random colors (reference random_colors)."""
    brightness = 1.0 if bright else 0.7
    hsv = [(i / N, 1, brightness) for i in range(N)]
    colors = list(map(lambda c: colorsys.hsv_to_rgb(*c), hsv))
    random.shuffle(colors)
    return colors


def apply_mask(image, mask, color, alpha=0.5):
    """Alpha-blend a binary mask onto an image (reference apply_mask)."""
    for c in range(3):
        image[:, :, c] = np.where(
            mask == 1,
            image[:, :, c] * (1 - alpha) + alpha * color[c] * 255,
            image[:, :, c])
    return image


def display_images(images, titles=None, cols=4, cmap=None, norm=None,
                   interpolation=None, save_path=None):
    plt = _plt()
    titles = titles if titles is not None else [""] * len(images)
    rows = len(images) // cols + 1
    plt.figure(figsize=(14, 14 * rows // cols))
    i = 1
    for image, title in zip(images, titles):
        plt.subplot(rows, cols, i)
        plt.title(title, fontsize=9)
        plt.axis('off')
        plt.imshow(image.astype(np.uint8), cmap=cmap, norm=norm,
                   interpolation=interpolation)
        i += 1
    if save_path:
        plt.savefig(save_path)
        plt.close()
    else:
        plt.show()


def display_instances(image, boxes, masks, class_ids, class_names,
                      scores=None, title="", figsize=(16, 16), ax=None,
                      show_mask=True, show_bbox=True, colors=None,
                      captions=None, save_path=None):
    """Draw boxes + masks + captions (reference display_instances).

    boxes: [N, (y1,x1,y2,x2)] pixel; masks: [H, W, N]; class_ids: [N].
    """
    plt = _plt()
    from matplotlib import patches

    N = boxes.shape[0]
    if not N:
        print("\n*** No instances to display *** \n")
    else:
        assert boxes.shape[0] == masks.shape[-1] == class_ids.shape[0]

    auto_show = False
    if not ax:
        _, ax = plt.subplots(1, figsize=figsize)
        auto_show = True

    colors = colors or random_colors(N)
    height, width = image.shape[:2]
    ax.set_ylim(height + 10, -10)
    ax.set_xlim(-10, width + 10)
    ax.axis('off')
    ax.set_title(title)

    masked_image = image.astype(np.uint32).copy()
    for i in range(N):
        color = colors[i]
        if not np.any(boxes[i]):
            continue
        y1, x1, y2, x2 = boxes[i]
        if show_bbox:
            p = patches.Rectangle((x1, y1), x2 - x1, y2 - y1, linewidth=2,
                                  alpha=0.7, linestyle="dashed",
                                  edgecolor=color, facecolor='none')
            ax.add_patch(p)

        if captions is None:
            class_id = class_ids[i]
            score = scores[i] if scores is not None else None
            label = class_names[class_id]
            caption = f"{label} {score:.3f}" if score else label
        else:
            caption = captions[i]
        ax.text(x1, y1 + 8, caption, color='w', size=11,
                backgroundcolor="none")

        mask = masks[:, :, i]
        if show_mask:
            masked_image = apply_mask(masked_image, mask, color)
        # Mask polygon outline.
        padded_mask = np.zeros(
            (mask.shape[0] + 2, mask.shape[1] + 2), dtype=np.uint8)
        padded_mask[1:-1, 1:-1] = mask
        for verts in find_contours(padded_mask, 0.5):
            verts = np.fliplr(verts) - 1
            p = patches.Polygon(verts, facecolor="none", edgecolor=color)
            ax.add_patch(p)
    ax.imshow(masked_image.astype(np.uint8))
    if save_path:
        plt.savefig(save_path, bbox_inches="tight")
        plt.close()
    elif auto_show:
        plt.show()
    return ax


def display_differences(image, gt_box, gt_class_id, gt_mask, pred_box,
                        pred_class_id, pred_score, pred_mask, class_names,
                        title="", ax=None, show_mask=True, show_box=True,
                        iou_threshold=0.5, score_threshold=0.5,
                        save_path=None):
    """GT (green) vs predictions (red/captioned) — reference parity."""
    gt_match, pred_match, overlaps = compute_matches(
        gt_box, gt_class_id, gt_mask, pred_box, pred_class_id, pred_score,
        pred_mask, iou_threshold=iou_threshold,
        score_threshold=score_threshold)
    colors = [(0, 1, 0, .8)] * len(gt_box) \
        + [(1, 0, 0, 1)] * len(pred_box)
    class_ids = np.concatenate([gt_class_id, pred_class_id])
    scores = np.concatenate([np.zeros([len(gt_box)]), pred_score])
    boxes = np.concatenate([gt_box, pred_box])
    masks = np.concatenate([gt_mask, pred_mask], axis=-1)
    captions = ["" for _ in range(len(gt_box))] + [
        "{:.2f} / {:.2f}".format(
            pred_score[i],
            (overlaps[i, int(pred_match[i])]
             if pred_match[i] > -1 else overlaps[i].max()))
        for i in range(len(pred_box))]
    title = title or "Ground Truth and Detections\n GT=green, pred=red, captions: score/IoU"
    return display_instances(
        image, boxes, masks, class_ids, class_names, scores, ax=ax,
        show_bbox=show_box, show_mask=show_mask, colors=colors,
        captions=captions, title=title, save_path=save_path)


def draw_rois(image, rois, refined_rois, mask, class_ids, class_names,
              limit=10, save_path=None):
    """Anchors/refined-roi debugging plot (reference draw_rois)."""
    plt = _plt()
    from matplotlib import patches

    masked_image = image.copy()
    ids = np.arange(rois.shape[0], dtype=np.int32)
    ids = np.random.choice(ids, limit,
                           replace=False) if ids.shape[0] > limit else ids
    fig, ax = plt.subplots(1, figsize=(12, 12))
    ax.set_ylim(image.shape[0] + 20, -20)
    ax.set_xlim(-50, image.shape[1] + 20)
    ax.axis('off')
    for i, id_ in enumerate(ids):
        color = np.random.rand(3)
        class_id = class_ids[id_]
        y1, x1, y2, x2 = rois[id_]
        p = patches.Rectangle((x1, y1), x2 - x1, y2 - y1, linewidth=2,
                              edgecolor=color if class_id else "gray",
                              facecolor='none', linestyle="dashed")
        ax.add_patch(p)
        if class_id:
            ry1, rx1, ry2, rx2 = refined_rois[id_]
            p = patches.Rectangle((rx1, ry1), rx2 - rx1, ry2 - ry1,
                                  linewidth=2, edgecolor=color,
                                  facecolor='none')
            ax.add_patch(p)
            ax.add_line(plt.matplotlib.lines.Line2D(
                [x1, rx1], [y1, ry1], color=color))
            label = class_names[class_id]
            ax.text(rx1, ry1 + 8, label, color='w', size=11,
                    backgroundcolor="none")
    ax.imshow(masked_image.astype(np.uint8))
    if save_path:
        plt.savefig(save_path)
        plt.close()


def draw_box(image, box, color):
    """In-place 2px box painting on a numpy image (reference draw_box)."""
    y1, x1, y2, x2 = box
    image[y1:y1 + 2, x1:x2] = color
    image[y2:y2 + 2, x1:x2] = color
    image[y1:y2, x1:x1 + 2] = color
    image[y1:y2, x2:x2 + 2] = color
    return image


def display_top_masks(image, mask, class_ids, class_names, limit=4,
                      save_path=None):
    """Largest masks per class (reference display_top_masks)."""
    to_display = []
    titles = []
    to_display.append(image)
    titles.append(f"H x W={image.shape[0]} x {image.shape[1]}")
    unique_class_ids = np.unique(class_ids)
    mask_area = [np.sum(mask[:, :, np.where(class_ids == i)[0]])
                 for i in unique_class_ids]
    top_ids = [v[0] for v in sorted(zip(unique_class_ids, mask_area),
                                    key=lambda r: r[1], reverse=True)
               if v[1] > 0]
    for i in range(limit):
        class_id = top_ids[i] if i < len(top_ids) else -1
        m = mask[:, :, np.where(class_ids == class_id)[0]]
        m = np.sum(m * np.arange(1, m.shape[-1] + 1), -1)
        to_display.append(m)
        titles.append(class_names[class_id] if class_id != -1 else "-")
    display_images(to_display, titles=titles, cols=limit + 1,
                   cmap="Blues_r", save_path=save_path)


def plot_precision_recall(AP, precisions, recalls, save_path=None):
    plt = _plt()
    _, ax = plt.subplots(1)
    ax.set_title(f"Precision-Recall Curve. AP@50 = {AP:.3f}")
    ax.set_ylim(0, 1.1)
    ax.set_xlim(0, 1.1)
    ax.plot(recalls, precisions)
    if save_path:
        plt.savefig(save_path)
        plt.close()


def plot_overlaps(gt_class_ids, pred_class_ids, pred_scores, overlaps,
                  class_names, threshold=0.5, save_path=None):
    """Grid of IoU overlaps pred x gt (reference plot_overlaps)."""
    plt = _plt()
    plt.figure(figsize=(12, 10))
    plt.imshow(overlaps, interpolation='nearest', cmap=plt.cm.Blues)
    plt.yticks(np.arange(len(pred_class_ids)),
               [f"{class_names[int(i)]} ({s:.2f})"
                for i, s in zip(pred_class_ids, pred_scores)])
    plt.xticks(np.arange(len(gt_class_ids)),
               [class_names[int(i)] for i in gt_class_ids], rotation=90)
    for i in range(overlaps.shape[0]):
        for j in range(overlaps.shape[1]):
            text = ""
            if overlaps[i, j] > threshold:
                text = "match" if gt_class_ids[j] == pred_class_ids[i] \
                    else "wrong"
            plt.text(j, i, f"{overlaps[i, j]:.3f}\n{text}",
                     horizontalalignment="center",
                     verticalalignment="center",
                     fontsize=9)
    plt.xlabel("Ground Truth")
    plt.ylabel("Predictions")
    if save_path:
        plt.savefig(save_path)
        plt.close()


def draw_boxes(image, boxes=None, refined_boxes=None, masks=None,
               captions=None, visibilities=None, title="", ax=None,
               save_path=None):
    """General-purpose box/mask drawing (reference draw_boxes)."""
    plt = _plt()
    from matplotlib import patches

    N = boxes.shape[0] if boxes is not None else \
        (refined_boxes.shape[0] if refined_boxes is not None else 0)
    auto_show = False
    if not ax:
        _, ax = plt.subplots(1, figsize=(12, 12))
        auto_show = True
    colors = random_colors(N)
    margin = image.shape[0] // 10
    ax.set_ylim(image.shape[0] + margin, -margin)
    ax.set_xlim(-margin, image.shape[1] + margin)
    ax.axis('off')
    ax.set_title(title)
    masked_image = image.astype(np.uint32).copy()
    for i in range(N):
        visibility = visibilities[i] if visibilities is not None else 1
        if visibility == 0:
            color, style, alpha = "gray", "dotted", 0.5
        elif visibility == 1:
            color, style, alpha = colors[i], "dotted", 1
        else:
            color, style, alpha = colors[i], "solid", 1
        if boxes is not None:
            if not np.any(boxes[i]):
                continue
            y1, x1, y2, x2 = boxes[i]
            p = patches.Rectangle((x1, y1), x2 - x1, y2 - y1, linewidth=2,
                                  alpha=alpha, linestyle=style,
                                  edgecolor=color, facecolor='none')
            ax.add_patch(p)
        if refined_boxes is not None and visibility > 0:
            ry1, rx1, ry2, rx2 = refined_boxes[i].astype(np.int32)
            p = patches.Rectangle((rx1, ry1), rx2 - rx1, ry2 - ry1,
                                  linewidth=2, edgecolor=color,
                                  facecolor='none')
            ax.add_patch(p)
            if boxes is not None:
                ax.add_line(plt.matplotlib.lines.Line2D(
                    [x1, rx1], [y1, ry1], color=color))
        if captions is not None:
            x = boxes[i][1] if boxes is not None else refined_boxes[i][1]
            y = boxes[i][0] if boxes is not None else refined_boxes[i][0]
            ax.text(x, y, captions[i], size=11, verticalalignment='top',
                    color='w', backgroundcolor="none")
        if masks is not None:
            mask = masks[:, :, i]
            masked_image = apply_mask(masked_image, mask, color)
    ax.imshow(masked_image.astype(np.uint8))
    if save_path:
        plt.savefig(save_path)
        plt.close()
    elif auto_show:
        plt.show()


def display_table(table):
    """Print a list-of-lists as a text table (reference display_table
    used IPython HTML; terminal-native here)."""
    widths = [max(len(str(row[i])) for row in table)
              for i in range(len(table[0]))]
    for row in table:
        print("  ".join(str(v).ljust(w) for v, w in zip(row, widths)))


def display_weight_stats(model):
    """Weight shape/min/max/std table (reference display_weight_stats).

    Accepts the MaskRCNN facade (reference signature) or a raw
    nn.Module."""
    net = getattr(model, "model", model)
    table = [["WEIGHT NAME", "SHAPE", "MIN", "MAX", "STD"]]
    for name, p in net.named_parameters():
        weight = p.detach().cpu().numpy()
        alert = ""
        if weight.min() == weight.max():
            alert = "*** dead?"
        if np.abs(weight).max() > 1000:
            alert += " *** overflow?"
        table.append([
            name + alert, str(tuple(weight.shape)),
            f"{weight.min():+9.4f}", f"{weight.max():+10.4f}",
            f"{weight.std():+9.4f}",
        ])
    display_table(table)
    return table
