"""URL resolution (parity: reference tests/test_fs_utils.py)."""
import pytest

from petastorm_amd.fs_utils import (get_filesystem_and_path_or_paths,
                                    normalize_dataset_url_or_urls,
                                    normalize_dir_url)


def test_normalize_dir_url():
    assert normalize_dir_url('file:///a/b/') == 'file:///a/b'
    with pytest.raises(ValueError):
        normalize_dir_url(123)


def test_normalize_url_list():
    assert normalize_dataset_url_or_urls(['file:///a/', 'file:///b']) == \
        ['file:///a', 'file:///b']
    with pytest.raises(ValueError):
        normalize_dataset_url_or_urls([])


def test_file_url_resolution(tmp_path):
    fs, path = get_filesystem_and_path_or_paths('file://' + str(tmp_path))
    assert path == str(tmp_path)
    assert fs.exists(str(tmp_path))


def test_bare_path_resolution(tmp_path):
    fs, path = get_filesystem_and_path_or_paths(str(tmp_path))
    assert path == str(tmp_path)


def test_mixed_schemes_raise():
    with pytest.raises(ValueError):
        get_filesystem_and_path_or_paths(['file:///a', 's3://bucket/b'])


def test_url_list_resolution(tmp_path):
    fs, paths = get_filesystem_and_path_or_paths(
        ['file://' + str(tmp_path), 'file://' + str(tmp_path)])
    assert paths == [str(tmp_path)] * 2
