"""Terminal plot of a logs.tsv column over steps.

Capability parity with the reference's examples/plot.py (gnuplot-style
terminal plots of training logs).

Usage: python examples/plot.py path/to/logs.tsv [column] [--width N] [--height N]
"""
import argparse
import csv


def load(path, column):
    xs, ys = [], []
    with open(path) as f:
        for row in csv.DictReader(f, delimiter="\t"):
            try:
                x = float(row.get("step", len(xs)))
                y = float(row[column])
            except (TypeError, ValueError, KeyError):
                continue
            xs.append(x)
            ys.append(y)
    return xs, ys


def plot(xs, ys, width=100, height=24, title=""):
    if not xs:
        print("(no data)")
        return
    xmin, xmax = min(xs), max(xs)
    ymin, ymax = min(ys), max(ys)
    if xmax == xmin:
        xmax += 1
    if ymax == ymin:
        ymax += 1
    grid = [[" "] * width for _ in range(height)]
    for x, y in zip(xs, ys):
        col = int((x - xmin) / (xmax - xmin) * (width - 1))
        row = int((y - ymin) / (ymax - ymin) * (height - 1))
        grid[height - 1 - row][col] = "*"
    print(title)
    for i, line in enumerate(grid):
        label = ymax - (ymax - ymin) * i / (height - 1)
        print("%12.2f |%s" % (label, "".join(line)))
    print(" " * 13 + "+" + "-" * width)
    print(" " * 14 + "%-.0f%s%.0f" % (xmin, " " * (width - 20), xmax))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("path")
    ap.add_argument("column", nargs="?", default="mean_episode_return")
    ap.add_argument("--width", type=int, default=100)
    ap.add_argument("--height", type=int, default=24)
    args = ap.parse_args()
    xs, ys = load(args.path, args.column)
    plot(xs, ys, args.width, args.height, title="%s (%s)" % (args.column, args.path))


if __name__ == "__main__":
    main()
