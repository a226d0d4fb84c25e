"""Convert any PIL-readable image to binary PPM (P6) for helmet_infer.

python tools/cpp_infer/to_ppm.py input.jpg [output.ppm]
"""

import sys

from PIL import Image

if __name__ == '__main__':
    src = sys.argv[1]
    dst = sys.argv[2] if len(sys.argv) > 2 else src.rsplit('.', 1)[0] + '.ppm'
    Image.open(src).convert('RGB').save(dst, format='PPM')
    print(dst)
