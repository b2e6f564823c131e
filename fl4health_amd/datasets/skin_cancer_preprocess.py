"""Raw-metadata preprocessor for the skin-cancer federation (capability of
reference fl4health/datasets/skin_cancer/preprocess_skin.py:1-332): turns each
public dataset's metadata CSV into a unified per-site JSON manifest whose rows
carry an image path plus one-hot labels in both the site's ORIGINAL diagnosis
space and the shared OFFICIAL 8-class space (MEL/NV/BCC/AK/BKL/DF/VASC/SCC).

Design differs from the reference (four near-identical per-dataset functions):
each site is a declarative ``SiteSpec`` — metadata location, image-path
template, diagnosis column, label map, native column list — and one engine
(`preprocess_site`) does the work. ISIC-2019 rows are already one-hot in the
CSV, so its spec uses `onehot_columns` instead of a label map; the HAM10000
CSV splits into two sites (rosendahl / vienna) on its `dataset` column.

Manifests feed `skin_cancer.load_skin_cancer_data` when real data is present.
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Callable

import pandas as pd

OFFICIAL_COLUMNS = ["MEL", "NV", "BCC", "AK", "BKL", "DF", "VASC", "SCC"]

HAM10000_LABEL_MAP = {
    "akiec": "AK",
    "bcc": "BCC",
    "bkl": "BKL",
    "df": "DF",
    "mel": "MEL",
    "nv": "NV",
    "vasc": "VASC",
}

PAD_UFES_20_LABEL_MAP = {
    "ACK": "AK",
    "BCC": "BCC",
    "MEL": "MEL",
    "NEV": "NV",
    "SCC": "SCC",
    "SEK": "BKL",
}

DERM7PT_LABEL_MAP = {
    "basal cell carcinoma": "BCC",
    "blue nevus": "NV",
    "clark nevus": "NV",
    "combined nevus": "NV",
    "congenital nevus": "NV",
    "dermal nevus": "NV",
    "dermatofibroma": "DF",
    "lentigo": "MISC",
    "melanoma": "MEL",
    "melanoma (0.76 to 1.5 mm)": "MEL",
    "melanoma (in situ)": "MEL",
    "melanoma (less than 0.76 mm)": "MEL",
    "melanoma (more than 1.5 mm)": "MEL",
    "melanoma metastasis": "MEL",
    "melanosis": "MISC",
    "miscellaneous": "MISC",
    "recurrent nevus": "NV",
    "reed or spitz nevus": "NV",
    "seborrheic keratosis": "BKL",
    "vascular lesion": "VASC",
}


@dataclass
class SiteSpec:
    """Everything the manifest engine needs to know about one federated site."""

    name: str  # manifest stem, e.g. "HAM_rosendahl"
    metadata: str  # CSV path relative to the dataset root
    image_dir: str  # joined with the image column to form img_path
    image_column: str  # CSV column holding the image id / filename
    image_suffix: str = ""  # appended to the image column value (".jpg" ...)
    diagnosis_column: str | None = None  # column mapped through label_map
    label_map: dict[str, str] = field(default_factory=dict)
    native_columns: list[str] = field(default_factory=list)
    onehot_columns: list[str] | None = None  # CSV already one-hot (ISIC-2019)
    row_filter: Callable[[pd.DataFrame], pd.DataFrame] | None = None


def save_to_json(data: dict[str, Any], path: str | Path) -> None:
    with open(path, "w", encoding="utf-8") as f:
        json.dump(data, f, indent="\t")


def one_hot(label: str, columns: list[str]) -> list[int]:
    """One-hot `label` against `columns`; unknown labels (e.g. MISC lesions
    outside the official 8) one-hot to all-zeros in that space."""
    return [1 if c == label else 0 for c in columns]


def preprocess_site(
    spec: SiteSpec, data_root: str | Path, official_columns: list[str] | None = None
) -> Path:
    """Build `<data_root>/<site>.json`: rows of {img_path, origin_labels
    (site-native one-hot), extended_labels (official-space one-hot)}."""
    official = official_columns or OFFICIAL_COLUMNS
    root = Path(data_root)
    df = pd.read_csv(root / spec.metadata)
    if spec.row_filter is not None:
        df = spec.row_filter(df)
    manifest: dict[str, Any] = {
        "columns": official,
        "original_columns": spec.onehot_columns or spec.native_columns,
        "data": [],
    }
    for _, row in df.iterrows():
        img = os.path.join(str(root), spec.image_dir, str(row[spec.image_column]) + spec.image_suffix)
        if spec.onehot_columns is not None:
            origin = [int(row[c]) for c in spec.onehot_columns]
            extended = [int(row[c]) if c in spec.onehot_columns else 0 for c in official]
        else:
            label = spec.label_map[row[spec.diagnosis_column]]
            origin = one_hot(label, spec.native_columns)
            extended = one_hot(label, official)
        manifest["data"].append(
            {"img_path": img, "origin_labels": origin, "extended_labels": extended}
        )
    out = root / f"{spec.name}.json"
    save_to_json(manifest, out)
    return out


def extract_isic_barcelona(data_root: str | Path) -> Path:
    """ISIC-2019's federated site is the Barcelona (BCN lesion-id) subset.
    Select it via the companion metadata CSV and write `ISIC_2019_core.csv`,
    which the ISIC SiteSpec then consumes as its metadata."""
    root = Path(data_root)
    gt = pd.read_csv(root / "ISIC_2019/ISIC_2019_Training_GroundTruth.csv")
    meta = pd.read_csv(root / "ISIC_2019/ISIC_2019_Training_Metadata.csv")
    bcn_ids = meta[meta["lesion_id"].fillna("").str.contains("BCN")]["image"]
    core = gt[gt["image"].isin(bcn_ids)].reset_index(drop=True)
    out = root / "ISIC_2019/ISIC_2019_core.csv"
    core.to_csv(out, index=False)
    return out


def default_site_specs() -> dict[str, SiteSpec]:
    """The four public datasets as six federated sites (ISIC counts once here;
    HAM10000 splits in two; reference preprocess_skin.py __main__:320-332)."""
    return {
        "ISIC_19_Barcelona": SiteSpec(
            name="ISIC_19_Barcelona",
            metadata="ISIC_2019/ISIC_2019_core.csv",
            image_dir="ISIC_2019/ISIC_2019_Training_Input",
            image_column="image",
            image_suffix=".jpg",
            onehot_columns=OFFICIAL_COLUMNS,
        ),
        "HAM_rosendahl": SiteSpec(
            name="HAM_rosendahl",
            metadata="HAM10000/HAM10000_metadata",
            image_dir="HAM10000",
            image_column="image_id",
            image_suffix=".jpg",
            diagnosis_column="dx",
            label_map=HAM10000_LABEL_MAP,
            native_columns=["MEL", "NV", "BCC", "AK", "BKL", "DF", "VASC"],
            row_filter=lambda df: df[df["dataset"] == "rosendahl"].reset_index(drop=True),
        ),
        "HAM_vienna": SiteSpec(
            name="HAM_vienna",
            metadata="HAM10000/HAM10000_metadata",
            image_dir="HAM10000",
            image_column="image_id",
            image_suffix=".jpg",
            diagnosis_column="dx",
            label_map=HAM10000_LABEL_MAP,
            native_columns=["MEL", "NV", "BCC", "AK", "BKL", "DF", "VASC"],
            row_filter=lambda df: df[df["dataset"] != "rosendahl"].reset_index(drop=True),
        ),
        "PAD_UFES_20": SiteSpec(
            name="PAD_UFES_20",
            metadata="PAD-UFES-20/metadata.csv",
            image_dir="PAD-UFES-20",
            image_column="img_id",
            diagnosis_column="diagnostic",
            label_map=PAD_UFES_20_LABEL_MAP,
            native_columns=["MEL", "NV", "BCC", "AK", "BKL", "SCC"],
        ),
        "Derm7pt": SiteSpec(
            name="Derm7pt",
            metadata="Derm7pt/meta/meta_core.csv",
            image_dir="Derm7pt/images",
            image_column="derm",
            diagnosis_column="diagnosis",
            label_map=DERM7PT_LABEL_MAP,
            native_columns=["MEL", "NV", "BCC", "BKL", "DF", "VASC"],
        ),
    }


def preprocess_all(data_root: str | Path, sites: list[str] | None = None) -> list[Path]:
    """Preprocess every requested site whose metadata exists under
    `data_root`; returns the manifests written. Missing sites are skipped
    (offline image: there is no downloaded data here)."""
    specs = default_site_specs()
    root = Path(data_root)
    written: list[Path] = []
    for name in sites or list(specs):
        spec = specs[name]
        if name == "ISIC_19_Barcelona" and (root / "ISIC_2019/ISIC_2019_Training_GroundTruth.csv").exists():
            extract_isic_barcelona(root)
        if not (root / spec.metadata).exists():
            continue
        written.append(preprocess_site(spec, root))
    return written
