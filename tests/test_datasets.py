

# ---------------------------------------------------------------------------
# skin-cancer raw-metadata preprocessor (reference preprocess_skin.py parity)
# ---------------------------------------------------------------------------

def _write_skin_fixture(root):
    """Synthetic metadata trees for all four datasets (six sites)."""
    import pandas as pd

    (root / "ISIC_2019").mkdir(parents=True)
    pd.DataFrame(
        {
            "image": ["ISIC_001", "ISIC_002", "ISIC_003"],
            "MEL": [1, 0, 0], "NV": [0, 1, 0], "BCC": [0, 0, 1],
            "AK": [0, 0, 0], "BKL": [0, 0, 0], "DF": [0, 0, 0],
            "VASC": [0, 0, 0], "SCC": [0, 0, 0], "UNK": [0, 0, 0],
        }
    ).to_csv(root / "ISIC_2019/ISIC_2019_Training_GroundTruth.csv", index=False)
    pd.DataFrame(
        {"image": ["ISIC_001", "ISIC_002", "ISIC_003"],
         "lesion_id": ["BCN_1", None, "HAM_9"]}
    ).to_csv(root / "ISIC_2019/ISIC_2019_Training_Metadata.csv", index=False)
    (root / "HAM10000").mkdir()
    pd.DataFrame(
        {"image_id": ["h1", "h2", "h3"], "dx": ["mel", "bkl", "akiec"],
         "dataset": ["rosendahl", "vienna", "vidir_modern"]}
    ).to_csv(root / "HAM10000/HAM10000_metadata", index=False)
    (root / "PAD-UFES-20").mkdir()
    pd.DataFrame(
        {"img_id": ["p1.png", "p2.png"], "diagnostic": ["ACK", "SEK"]}
    ).to_csv(root / "PAD-UFES-20/metadata.csv", index=False)
    (root / "Derm7pt/meta").mkdir(parents=True)
    pd.DataFrame(
        {"derm": ["d1.jpg", "d2.jpg", "d3.jpg"],
         "diagnosis": ["melanoma (in situ)", "clark nevus", "lentigo"]}
    ).to_csv(root / "Derm7pt/meta/meta_core.csv", index=False)


def test_skin_cancer_preprocess_all_sites(tmp_path):
    import json

    from fl4health_amd.datasets.skin_cancer_preprocess import (
        OFFICIAL_COLUMNS,
        preprocess_all,
    )

    _write_skin_fixture(tmp_path)
    manifests = {p.stem: p for p in preprocess_all(tmp_path)}
    assert set(manifests) == {
        "ISIC_19_Barcelona", "HAM_rosendahl", "HAM_vienna", "PAD_UFES_20", "Derm7pt"
    }
    # ISIC: only the BCN lesion survives the Barcelona filter, one-hot intact
    isic = json.loads(manifests["ISIC_19_Barcelona"].read_text())
    assert len(isic["data"]) == 1
    assert isic["data"][0]["img_path"].endswith("ISIC_001.jpg")
    assert isic["data"][0]["extended_labels"] == [1, 0, 0, 0, 0, 0, 0, 0]
    # HAM splits on the dataset column: rosendahl=1 row, vienna=2 rows
    ros = json.loads(manifests["HAM_rosendahl"].read_text())
    vie = json.loads(manifests["HAM_vienna"].read_text())
    assert len(ros["data"]) == 1 and len(vie["data"]) == 2
    assert ros["data"][0]["extended_labels"][OFFICIAL_COLUMNS.index("MEL")] == 1
    # label remap: HAM akiec->AK; native space is 7-wide, official 8-wide
    akiec = vie["data"][1]
    assert akiec["extended_labels"][OFFICIAL_COLUMNS.index("AK")] == 1
    assert len(akiec["origin_labels"]) == 7 and len(akiec["extended_labels"]) == 8
    # PAD: ACK->AK, SEK->BKL
    pad = json.loads(manifests["PAD_UFES_20"].read_text())
    assert pad["data"][0]["extended_labels"][OFFICIAL_COLUMNS.index("AK")] == 1
    assert pad["data"][1]["extended_labels"][OFFICIAL_COLUMNS.index("BKL")] == 1
    assert pad["data"][0]["img_path"].endswith("p1.png")  # no suffix appended
    # Derm7pt: melanoma variants->MEL, nevus variants->NV, MISC -> all-zero
    derm = json.loads(manifests["Derm7pt"].read_text())
    assert derm["data"][0]["extended_labels"][OFFICIAL_COLUMNS.index("MEL")] == 1
    assert derm["data"][1]["extended_labels"][OFFICIAL_COLUMNS.index("NV")] == 1
    assert sum(derm["data"][2]["extended_labels"]) == 0  # lentigo = MISC


def test_skin_cancer_preprocess_skips_missing_sites(tmp_path):
    from fl4health_amd.datasets.skin_cancer_preprocess import preprocess_all

    assert preprocess_all(tmp_path) == []  # empty root: nothing written, no raise
