"""Amino-acid constants for the all-atom geometry.

Standard public biochemistry data (IUPAC atom nomenclature, side-chain
chi-angle definitions) in the AlphaFold atom37 convention — covers the
tables the torsion/frames code needs from the reference's
residue_constants.py (ppfleetx/models/protein_folding/
residue_constants.py, itself standard AlphaFold data).
"""

from __future__ import annotations

# 20 restypes, alphabetical by 3-letter code (the AlphaFold order)
restypes = [
    "A", "R", "N", "D", "C", "Q", "E", "G", "H", "I",
    "L", "K", "M", "F", "P", "S", "T", "W", "Y", "V",
]
restype_order = {r: i for i, r in enumerate(restypes)}
restype_num = len(restypes)  # 20; index 20 = UNK

restype_1to3 = {
    "A": "ALA", "R": "ARG", "N": "ASN", "D": "ASP", "C": "CYS",
    "Q": "GLN", "E": "GLU", "G": "GLY", "H": "HIS", "I": "ILE",
    "L": "LEU", "K": "LYS", "M": "MET", "F": "PHE", "P": "PRO",
    "S": "SER", "T": "THR", "W": "TRP", "Y": "TYR", "V": "VAL",
}
restype_3to1 = {v: k for k, v in restype_1to3.items()}

# the 37 heavy-atom types (atom37 convention), fixed order
atom_types = [
    "N", "CA", "C", "CB", "O", "CG", "CG1", "CG2", "OG", "OG1", "SG",
    "CD", "CD1", "CD2", "ND1", "ND2", "OD1", "OD2", "SD", "CE", "CE1",
    "CE2", "CE3", "NE", "NE1", "NE2", "OE1", "OE2", "CH2", "NH1", "NH2",
    "OH", "CZ", "CZ2", "CZ3", "NZ", "OXT",
]
atom_order = {a: i for i, a in enumerate(atom_types)}
atom_type_num = len(atom_types)  # 37

# side-chain chi angles: 4 defining atoms each, per 3-letter restype
chi_angles_atoms = {
    "ALA": [],
    "ARG": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD"],
            ["CB", "CG", "CD", "NE"], ["CG", "CD", "NE", "CZ"]],
    "ASN": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "OD1"]],
    "ASP": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "OD1"]],
    "CYS": [["N", "CA", "CB", "SG"]],
    "GLN": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD"],
            ["CB", "CG", "CD", "OE1"]],
    "GLU": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD"],
            ["CB", "CG", "CD", "OE1"]],
    "GLY": [],
    "HIS": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "ND1"]],
    "ILE": [["N", "CA", "CB", "CG1"], ["CA", "CB", "CG1", "CD1"]],
    "LEU": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD1"]],
    "LYS": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD"],
            ["CB", "CG", "CD", "CE"], ["CG", "CD", "CE", "NZ"]],
    "MET": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "SD"],
            ["CB", "CG", "SD", "CE"]],
    "PHE": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD1"]],
    "PRO": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD"]],
    "SER": [["N", "CA", "CB", "OG"]],
    "THR": [["N", "CA", "CB", "OG1"]],
    "TRP": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD1"]],
    "TYR": [["N", "CA", "CB", "CG"], ["CA", "CB", "CG", "CD1"]],
    "VAL": [["N", "CA", "CB", "CG1"]],
}

# chi_angles_mask[restype][chi] = chi exists for that restype
chi_angles_mask = [
    [1.0 if c < len(chi_angles_atoms[restype_1to3[r]]) else 0.0
     for c in range(4)]
    for r in restypes
] + [[0.0, 0.0, 0.0, 0.0]]  # UNK

# chis that are pi-periodic (terminal group 180-degree symmetric):
# ASP chi2, GLU chi3, PHE chi2, TYR chi2
chi_pi_periodic = []
for r in restypes:
    r3name = restype_1to3[r]
    row = [0.0, 0.0, 0.0, 0.0]
    if r3name == "ASP":
        row[1] = 1.0
    elif r3name == "GLU":
        row[2] = 1.0
    elif r3name in ("PHE", "TYR"):
        row[1] = 1.0
    chi_pi_periodic.append(row)
chi_pi_periodic.append([0.0, 0.0, 0.0, 0.0])  # UNK

# per-restype atom37 existence mask
restype_name_to_atoms = {
    "ALA": ["N", "CA", "C", "CB", "O"],
    "ARG": ["N", "CA", "C", "CB", "O", "CG", "CD", "NE", "NH1", "NH2", "CZ"],
    "ASN": ["N", "CA", "C", "CB", "O", "CG", "ND2", "OD1"],
    "ASP": ["N", "CA", "C", "CB", "O", "CG", "OD1", "OD2"],
    "CYS": ["N", "CA", "C", "CB", "O", "SG"],
    "GLN": ["N", "CA", "C", "CB", "O", "CG", "CD", "NE2", "OE1"],
    "GLU": ["N", "CA", "C", "CB", "O", "CG", "CD", "OE1", "OE2"],
    "GLY": ["N", "CA", "C", "O"],
    "HIS": ["N", "CA", "C", "CB", "O", "CG", "CD2", "ND1", "CE1", "NE2"],
    "ILE": ["N", "CA", "C", "CB", "O", "CG1", "CG2", "CD1"],
    "LEU": ["N", "CA", "C", "CB", "O", "CG", "CD1", "CD2"],
    "LYS": ["N", "CA", "C", "CB", "O", "CG", "CD", "CE", "NZ"],
    "MET": ["N", "CA", "C", "CB", "O", "CG", "SD", "CE"],
    "PHE": ["N", "CA", "C", "CB", "O", "CG", "CD1", "CD2", "CE1", "CE2",
            "CZ"],
    "PRO": ["N", "CA", "C", "CB", "O", "CG", "CD"],
    "SER": ["N", "CA", "C", "CB", "O", "OG"],
    "THR": ["N", "CA", "C", "CB", "O", "CG2", "OG1"],
    "TRP": ["N", "CA", "C", "CB", "O", "CG", "CD1", "CD2", "CE2", "CE3",
            "NE1", "CH2", "CZ2", "CZ3"],
    "TYR": ["N", "CA", "C", "CB", "O", "CG", "CD1", "CD2", "CE1", "CE2",
            "OH", "CZ"],
    "VAL": ["N", "CA", "C", "CB", "O", "CG1", "CG2"],
}

restype_atom37_mask = [
    [1.0 if a in restype_name_to_atoms[restype_1to3[r]] else 0.0
     for a in atom_types]
    for r in restypes
] + [[0.0] * atom_type_num]  # UNK
