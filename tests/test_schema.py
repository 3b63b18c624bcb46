import pyarrow as pa
import pytest

from fugue_amd.schema import Schema, SchemaError, expression_to_schema, to_pa_datatype


def test_parse_simple():
    s = Schema("a:int,b:str")
    assert s.names == ["a", "b"]
    assert s.types == [pa.int32(), pa.string()]
    assert str(s) == "a:int,b:str"


def test_parse_all_types():
    s = Schema(
        "a:bool,b:byte,c:short,d:int,e:long,f:float,g:double,h:str,"
        "i:bytes,j:date,k:datetime,l:decimal(5,2)"
    )
    assert s["e"].type == pa.int64()
    assert s["k"].type == pa.timestamp("us")
    assert s["l"].type == pa.decimal128(5, 2)


def test_parse_nested():
    s = Schema("a:[int],b:{x:long,y:[str]},c:<str,int>")
    assert s["a"].type == pa.list_(pa.int32())
    assert s["b"].type == pa.struct([pa.field("x", pa.int64()), pa.field("y", pa.list_(pa.string()))])
    assert s["c"].type == pa.map_(pa.string(), pa.int32())
    # round trip
    assert Schema(str(s)) == s


def test_construct_variants():
    assert Schema(a=int, b=str) == "a:long,b:str"
    assert Schema([("a", "int"), ("b", pa.string())]) == "a:int,b:str"
    assert Schema(dict(a="int", b="str")) == "a:int,b:str"
    assert Schema("a:int", "b:str") == "a:int,b:str"
    assert Schema(Schema("a:int"), "b:str") == "a:int,b:str"


def test_errors():
    with pytest.raises(Exception):
        Schema("a:int,a:str")
    with pytest.raises(Exception):
        Schema("a:unknowntype")
    with pytest.raises(Exception):
        Schema(":int")


def test_ops():
    s = Schema("a:int,b:str,c:double")
    assert "a" in s
    assert "a:int" in s
    assert "a:str" not in s
    assert ["a", "b"] in s
    assert s.extract(["c", "a"]) == "c:double,a:int"
    assert s.exclude("b") == "a:int,c:double"
    assert s - "c:double" == "a:int,b:str"
    assert s + "d:bool" == "a:int,b:str,c:double,d:bool"
    assert s.rename({"a": "aa"}) == "aa:int,b:str,c:double"
    assert s.alter("a:long") == "a:long,b:str,c:double"
    assert s.union("b:str,d:int") == "a:int,b:str,c:double,d:int"
    assert s.intersect("c:double,b:str") == "b:str,c:double"


def test_uuid_stable():
    assert Schema("a:int").__uuid__() == Schema("a:int").__uuid__()
    assert Schema("a:int").__uuid__() != Schema("a:long").__uuid__()


def test_to_pa_datatype():
    assert to_pa_datatype("int") == pa.int32()
    assert to_pa_datatype(int) == pa.int64()
    assert to_pa_datatype("[long]") == pa.list_(pa.int64())
    assert to_pa_datatype(pa.int8()) == pa.int8()
