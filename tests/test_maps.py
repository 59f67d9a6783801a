"""Map type and functions (MapColumn offsets+keys+values layout)."""
import pytest

import sail_amd


@pytest.fixture()
def s():
    return sail_amd.SessionContext(device="cpu")


def test_map_construction(s):
    assert s.sql("SELECT map('a', 1, 'b', 2)").collect() == [({"a": 1, "b": 2},)]
    assert s.sql("SELECT map_from_arrays(array('x','y'), array(1,2))").collect() == [
        ({"x": 1, "y": 2},)]


def test_map_keys_values_size(s):
    rows = s.sql("SELECT map_keys(map('a',1)), map_values(map('a',1)), "
                 "size(map('a',1,'b',2))").collect()
    assert rows == [(["a"], [1], 2)]


def test_element_at_and_subscript(s):
    rows = s.sql("SELECT element_at(map('a',1,'b',2), 'b'), "
                 "element_at(map('a',1), 'z'), map('a',1)['a']").collect()
    assert rows == [(2, None, 1)]


def test_map_contains_key(s):
    rows = s.sql("SELECT map_contains_key(map('a',1), 'a'), "
                 "map_contains_key(map('a',1), 'z')").collect()
    assert rows == [(True, False)]


def test_per_row_key_lookup(s):
    s.create_dataframe({"k": ["p", "q"], "v": [1, 2]}, name="t")
    rows = s.sql("SELECT map(k, v)[k], map('p', 10, 'q', 20)[k] FROM t ORDER BY v").collect()
    assert rows == [(1, 10), (2, 20)]


def test_array_subscript_zero_based(s):
    assert s.sql("SELECT array(10, 20, 30)[1], array(10)[5]").collect() == [(20, None)]


def test_struct_roundtrip(s):
    s.create_dataframe({"a": [1, 2], "b": ["x", "y"]}, name="st")
    rows = s.sql("SELECT struct(a, b) FROM st ORDER BY a").collect()
    assert rows == [({"a": 1, "b": "x"},), ({"a": 2, "b": "y"},)]
    rows = s.sql("SELECT named_struct('p', a, 'q', b) FROM st ORDER BY a").collect()
    assert rows == [({"p": 1, "q": "x"},), ({"p": 2, "q": "y"},)]


def test_struct_field_access(s):
    s.create_dataframe({"a": [1, 2], "b": ["x", "y"]}, name="st2")
    assert s.sql("SELECT struct(a, b).b FROM st2 ORDER BY a").collect() == [("x",), ("y",)]
    assert s.sql("SELECT get_field(struct(a, b), 'a') + 10 FROM st2 ORDER BY a").collect() == [
        (11,), (12,)]
