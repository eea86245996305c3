#!/usr/bin/env python3
"""Annotation converters: voc<->yolo<->coco (reference: others/label_convert/ 8 scripts, folded into one CLI)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse
import json
import xml.etree.ElementTree as ET


def voc_to_yolo(xml_file, class_names):
    """VOC xml -> yolo txt lines (cls cx cy w h, normalized)."""
    tree = ET.parse(xml_file)
    size = tree.find("size")
    W = float(size.find("width").text)
    H = float(size.find("height").text)
    lines = []
    for obj in tree.findall("object"):
        name = obj.find("name").text
        if name not in class_names:
            continue
        cls = class_names.index(name)
        bb = obj.find("bndbox")
        x1, y1, x2, y2 = (float(bb.find(k).text)
                          for k in ("xmin", "ymin", "xmax", "ymax"))
        lines.append(f"{cls} {(x1+x2)/2/W:.6f} {(y1+y2)/2/H:.6f} "
                     f"{(x2-x1)/W:.6f} {(y2-y1)/H:.6f}")
    return lines


def yolo_to_voc_boxes(txt_lines, img_w, img_h):
    """yolo txt lines -> list of (cls, x1, y1, x2, y2) pixels."""
    out = []
    for line in txt_lines:
        c, cx, cy, w, h = (float(v) for v in line.split())
        out.append((int(c), (cx - w / 2) * img_w, (cy - h / 2) * img_h,
                    (cx + w / 2) * img_w, (cy + h / 2) * img_h))
    return out


def voc_to_coco(xml_files, class_names, out_json):
    """VOC xml set -> one COCO instances json."""
    images, annotations = [], []
    ann_id = 1
    for img_id, xf in enumerate(xml_files, 1):
        tree = ET.parse(xf)
        size = tree.find("size")
        W = int(size.find("width").text)
        H = int(size.find("height").text)
        fname = tree.find("filename").text
        images.append({"id": img_id, "file_name": fname,
                       "width": W, "height": H})
        for obj in tree.findall("object"):
            name = obj.find("name").text
            if name not in class_names:
                continue
            bb = obj.find("bndbox")
            x1, y1, x2, y2 = (float(bb.find(k).text)
                              for k in ("xmin", "ymin", "xmax", "ymax"))
            annotations.append({
                "id": ann_id, "image_id": img_id,
                "category_id": class_names.index(name) + 1,
                "bbox": [x1, y1, x2 - x1, y2 - y1],
                "area": (x2 - x1) * (y2 - y1), "iscrowd": 0})
            ann_id += 1
    coco = {"images": images, "annotations": annotations,
            "categories": [{"id": i + 1, "name": n}
                           for i, n in enumerate(class_names)]}
    with open(out_json, "w") as f:
        json.dump(coco, f)
    return coco


def yolo_to_voc_xml(txt_file, class_names, img_w, img_h, filename="img.jpg"):
    """yolo txt -> VOC annotation XML string."""
    from xml.dom import minidom
    from xml.etree.ElementTree import Element, SubElement, tostring

    root = Element("annotation")
    SubElement(root, "filename").text = filename
    size = SubElement(root, "size")
    SubElement(size, "width").text = str(img_w)
    SubElement(size, "height").text = str(img_h)
    SubElement(size, "depth").text = "3"
    with open(txt_file) as f:
        lines = [l for l in f if l.strip()]
    for cls, x1, y1, x2, y2 in yolo_to_voc_boxes(lines, img_w, img_h):
        obj = SubElement(root, "object")
        SubElement(obj, "name").text = class_names[cls]
        SubElement(obj, "difficult").text = "0"
        bb = SubElement(obj, "bndbox")
        SubElement(bb, "xmin").text = str(int(round(x1)))
        SubElement(bb, "ymin").text = str(int(round(y1)))
        SubElement(bb, "xmax").text = str(int(round(x2)))
        SubElement(bb, "ymax").text = str(int(round(y2)))
    return minidom.parseString(tostring(root)).toprettyxml(indent="  ")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("mode", choices=["voc2yolo", "voc2coco", "yolo2voc"])
    p.add_argument("inputs", nargs="+")
    p.add_argument("--classes", nargs="+", required=True)
    p.add_argument("--out", default="out.json")
    p.add_argument("--img-size", type=int, nargs=2, default=[640, 640],
                   help="width height (yolo2voc)")
    args = p.parse_args()
    if args.mode == "voc2yolo":
        for xf in args.inputs:
            print("\n".join(voc_to_yolo(xf, args.classes)))
    elif args.mode == "yolo2voc":
        for tf in args.inputs:
            print(yolo_to_voc_xml(tf, args.classes, *args.img_size))
    else:
        voc_to_coco(args.inputs, args.classes, args.out)
        print(f"wrote {args.out}")
